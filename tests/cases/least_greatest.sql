CREATE TABLE lg (ts TIMESTAMP TIME INDEX, host STRING PRIMARY KEY, a DOUBLE, b DOUBLE);
INSERT INTO lg VALUES (1000,'x',1,9),(2000,'x',7,3),(3000,'x',5,5);
SELECT ts, least(a,b), greatest(a,b) FROM lg ORDER BY ts;
SELECT greatest(least(a,b), 4) FROM lg ORDER BY ts;
