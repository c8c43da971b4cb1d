CREATE TABLE tzq (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO tzq (h, ts, v) VALUES ('a', 1451678400000, 1.0), ('a', 1451707200000, 2.0);
SELECT date_trunc('day', ts) AS d, count(*) AS c FROM tzq GROUP BY d ORDER BY d;
SET time_zone = '+08:00';
SELECT date_trunc('day', ts) AS d, count(*) AS c FROM tzq GROUP BY d ORDER BY d;
SHOW VARIABLES LIKE 'time_zone'
