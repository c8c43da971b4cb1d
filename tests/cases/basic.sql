CREATE TABLE g (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO g (h, ts, v) VALUES ('a', 1000, 1.0), ('b', 2000, 2.5), ('a', 3000, 3.0);
SELECT h, ts, v FROM g ORDER BY ts;
SELECT count(*), sum(v), min(v), max(v), avg(v) FROM g;
SELECT h, count(*) FROM g GROUP BY h ORDER BY h;
SELECT h, v FROM g WHERE v > 1.5 ORDER BY v;
SELECT h FROM g WHERE ts BETWEEN 1500 AND 2500;
