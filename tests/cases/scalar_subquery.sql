CREATE TABLE ssq (ts TIMESTAMP TIME INDEX, host STRING PRIMARY KEY, v DOUBLE);
INSERT INTO ssq VALUES (1000,'a',1),(2000,'b',5),(3000,'c',9);
SELECT host, v FROM ssq WHERE v > (SELECT avg(v) FROM ssq) ORDER BY host;
SELECT host, v - (SELECT min(v) FROM ssq) FROM ssq ORDER BY host;
