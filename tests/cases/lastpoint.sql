CREATE TABLE lp (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO lp (h, ts, v) VALUES ('a',1,1.0),('a',5,5.0),('a',3,3.0),('b',2,20.0),('b',9,90.0);
SELECT h, last_value(v) FROM lp GROUP BY h ORDER BY h;
SELECT h, max(ts) AS latest FROM lp GROUP BY h ORDER BY h
