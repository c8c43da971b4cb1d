CREATE TABLE er (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
SELECT count(*) FROM er;
SELECT h, v FROM er WHERE v > 100;
SELECT h, max(v) FROM er GROUP BY h;
INSERT INTO er (h, ts, v) VALUES ('a', 1, 1.0);
SELECT count(*) FROM er WHERE h = 'nope';
SELECT sum(v) FROM er WHERE h = 'nope'
