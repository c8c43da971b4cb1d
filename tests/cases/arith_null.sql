CREATE TABLE an (ts TIMESTAMP TIME INDEX, host STRING PRIMARY KEY, a DOUBLE, b DOUBLE);
INSERT INTO an VALUES (1000,'x',10,2),(2000,'x',5,0),(3000,'x',NULL,4);
SELECT ts, a + b, a - b, a * b FROM an ORDER BY ts;
SELECT ts, a / b FROM an ORDER BY ts;
SELECT sum(a), avg(b), count(a) FROM an;
