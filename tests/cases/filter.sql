CREATE TABLE f1 (h STRING, r STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h, r));
INSERT INTO f1 (h, r, ts, v) VALUES ('a','east',1,1.0),('b','west',2,2.0),('c','east',3,3.0),('d','west',4,4.0);
SELECT h FROM f1 WHERE v > 1.5 AND v < 3.5 ORDER BY h;
SELECT h FROM f1 WHERE r = 'east' ORDER BY h;
SELECT h FROM f1 WHERE r != 'east' ORDER BY h;
SELECT h FROM f1 WHERE h IN ('a', 'd') ORDER BY h;
SELECT h FROM f1 WHERE h NOT IN ('a', 'd') ORDER BY h;
SELECT h FROM f1 WHERE v BETWEEN 2 AND 3 ORDER BY h;
SELECT h FROM f1 WHERE h LIKE 'a%' OR h LIKE '%d' ORDER BY h;
SELECT h FROM f1 WHERE ts >= 2 AND ts <= 3 ORDER BY h
