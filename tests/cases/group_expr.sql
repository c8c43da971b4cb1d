CREATE TABLE ge (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO ge (h, ts, v) VALUES ('a',1000,1.0),('a',61000,2.0),('b',1000,3.0),('b',121000,4.0);
SELECT date_trunc('minute', ts) AS m, sum(v) AS s FROM ge GROUP BY m ORDER BY m;
SELECT h, date_trunc('minute', ts) AS m, max(v) AS mx FROM ge GROUP BY h, m ORDER BY h, m
