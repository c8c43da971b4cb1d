CREATE TABLE mp (ts TIMESTAMP TIME INDEX, dc STRING, host STRING, v DOUBLE, PRIMARY KEY (dc, host));
INSERT INTO mp VALUES (1000,'us','a',1),(2000,'us','b',2),(3000,'eu','a',3),(4000,'eu','c',4);
SELECT dc, host, v FROM mp ORDER BY dc, host;
SELECT dc, sum(v) FROM mp GROUP BY dc ORDER BY dc;
SELECT dc, host, max(v) FROM mp GROUP BY dc, host ORDER BY dc, host;
SELECT host, count(*) FROM mp WHERE dc = 'us' GROUP BY host ORDER BY host;
