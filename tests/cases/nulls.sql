CREATE TABLE n1 (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, w DOUBLE, PRIMARY KEY (h));
INSERT INTO n1 (h, ts, v, w) VALUES ('a',1,1.0,NULL),('b',2,NULL,2.0),('c',3,3.0,3.0);
SELECT h FROM n1 WHERE v IS NULL;
SELECT h FROM n1 WHERE v IS NOT NULL ORDER BY h;
SELECT count(v) AS cv, count(w) AS cw, count(*) AS call FROM n1;
SELECT sum(v) AS sv, avg(w) AS aw FROM n1;
SELECT h, coalesce(v, w, 0.0) AS c FROM n1 ORDER BY h
