CREATE TABLE hv (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h)) WITH ('append_mode'='true');
INSERT INTO hv (h, ts, v) VALUES ('a',1,1.0),('a',2,2.0),('b',3,10.0),('c',4,5.0);
SELECT h, sum(v) AS s FROM hv GROUP BY h HAVING sum(v) >= 5 ORDER BY h;
SELECT h, count(*) AS c FROM hv GROUP BY h HAVING count(*) > 1;
SELECT h, avg(v) AS a FROM hv GROUP BY h HAVING avg(v) < 2 AND count(*) = 2
