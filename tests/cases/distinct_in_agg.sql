CREATE TABLE da (h STRING, r STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h, r)) WITH ('append_mode'='true');
INSERT INTO da (h, r, ts, v) VALUES ('a','e',1,1.0),('a','w',2,1.0),('b','e',3,2.0),('b','e',4,3.0);
SELECT count(DISTINCT h) FROM da;
SELECT count(DISTINCT r) FROM da;
SELECT h, count(DISTINCT r) FROM da GROUP BY h ORDER BY h
