CREATE TABLE flv (ts TIMESTAMP TIME INDEX, host STRING PRIMARY KEY, v DOUBLE);
INSERT INTO flv VALUES (1000,'a',1),(3000,'a',3),(2000,'a',2),(1000,'b',10),(4000,'b',40);
SELECT host, last_value(v ORDER BY ts) FROM flv GROUP BY host ORDER BY host;
SELECT host, first_value(v ORDER BY ts) FROM flv GROUP BY host ORDER BY host;
