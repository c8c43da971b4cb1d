CREATE TABLE pq (ts TIMESTAMP TIME INDEX, dc STRING, h STRING, v DOUBLE, PRIMARY KEY (dc, h)) PARTITION ON COLUMNS (dc) (dc < 'g', dc >= 'g' AND dc < 'p', dc >= 'p');
INSERT INTO pq VALUES (1000,'ams','a',1),(2000,'lhr','b',2),(3000,'sfo','c',3),(4000,'ams','d',4);
SELECT dc, h, v FROM pq ORDER BY dc, h;
SELECT dc, sum(v) FROM pq GROUP BY dc ORDER BY dc;
SELECT count(*) FROM pq WHERE dc = 'ams';
SHOW CREATE TABLE pq;
