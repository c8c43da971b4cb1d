CREATE TABLE df (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO df VALUES (1000,'a',1),(61000,'a',2),(3601000,'a',3),(86401000,'a',4);
SELECT date_trunc('minute', ts), sum(v) FROM df GROUP BY date_trunc('minute', ts) ORDER BY 1;
SELECT date_trunc('hour', ts), count(*) FROM df GROUP BY date_trunc('hour', ts) ORDER BY 1;
SELECT date_bin(INTERVAL '1 hour', ts), max(v) FROM df GROUP BY date_bin(INTERVAL '1 hour', ts) ORDER BY 1;
