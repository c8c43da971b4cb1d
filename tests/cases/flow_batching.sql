CREATE TABLE fsrc (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
CREATE FLOW fgold SINK TO fsink AS SELECT h, date_bin('1 minute', ts) AS m, sum(v) AS s FROM fsrc GROUP BY h, m;
INSERT INTO fsrc (h, ts, v) VALUES ('a', 1000, 1.0), ('a', 2000, 2.0), ('b', 61000, 5.0);
ADMIN flow_tick();
SELECT h, m, s FROM fsink ORDER BY h
