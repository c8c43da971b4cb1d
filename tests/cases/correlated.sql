CREATE TABLE co (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h)) WITH ('append_mode'='true');
INSERT INTO co (h, ts, v) VALUES ('a',1,1.0),('a',2,3.0),('b',1,10.0),('b',2,20.0);
SELECT h, v FROM co t WHERE v > (SELECT avg(v) FROM co t2 WHERE t2.h = t.h) ORDER BY h;
SELECT count(*) FROM co t WHERE v = (SELECT max(v) FROM co t2 WHERE t2.h = t.h)
