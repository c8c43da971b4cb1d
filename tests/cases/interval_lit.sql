CREATE TABLE iv (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO iv (h, ts, v) VALUES ('a',0,1.0),('a',30000,2.0),('a',90000,3.0);
SELECT date_bin('1 minute', ts) AS m, count(*) AS c FROM iv GROUP BY m ORDER BY m;
SELECT date_bin('30 seconds', ts) AS m, sum(v) AS s FROM iv GROUP BY m ORDER BY m
