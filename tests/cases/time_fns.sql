CREATE TABLE tf (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO tf (h, ts, v) VALUES ('a', 1451606400000, 1.0), ('a', 1451610000000, 2.0), ('a', 1451692800000, 3.0);
SELECT date_trunc('hour', ts) AS hr, sum(v) AS s FROM tf GROUP BY hr ORDER BY hr;
SELECT date_trunc('day', ts) AS d, count(*) AS c FROM tf GROUP BY d ORDER BY d;
SELECT h, max(v) FROM tf WHERE ts >= 1451606400000 AND ts < 1451620000000 GROUP BY h
