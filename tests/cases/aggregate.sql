CREATE TABLE a1 (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO a1 (h, ts, v) VALUES ('x',1,1.0),('x',2,2.0),('y',1,10.0),('y',2,20.0),('y',3,NULL);
SELECT count(*) FROM a1;
SELECT count(v) FROM a1;
SELECT h, count(*) AS c, sum(v) AS s, avg(v) AS a, min(v) AS lo, max(v) AS hi FROM a1 GROUP BY h ORDER BY h;
SELECT sum(v) + count(*) AS mixed FROM a1;
SELECT h FROM a1 GROUP BY h HAVING sum(v) > 5 ORDER BY h;
SELECT count(DISTINCT h) FROM a1
