CREATE TABLE rw (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO rw (h, ts, v) VALUES ('a', 0, 1.0), ('a', 5000, 2.0), ('a', 10000, 4.0), ('a', 15000, 8.0), ('b', 0, 10.0), ('b', 10000, 20.0);
SELECT ts, h, sum(v) RANGE '10s' AS s FROM rw ALIGN '10s' ORDER BY h, ts;
SELECT ts, h, max(v) RANGE '20s' AS m FROM rw ALIGN '10s' ORDER BY h, ts;
SELECT ts, avg(v) RANGE '10s' AS a FROM rw ALIGN '10s' BY () ORDER BY ts;
SELECT ts, h, min(v) RANGE '5s' FILL PREV AS mn FROM rw WHERE h = 'b' ALIGN '5s' ORDER BY ts;
SELECT h, ts, row_number() OVER (PARTITION BY h ORDER BY ts) AS rn FROM rw ORDER BY h, ts;
SELECT h, ts, v - lag(v) OVER (PARTITION BY h ORDER BY ts) AS d FROM rw ORDER BY h, ts;
SELECT h, v / sum(v) OVER (PARTITION BY h) AS frac FROM rw ORDER BY h, v;
