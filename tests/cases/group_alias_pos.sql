CREATE TABLE gap (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO gap VALUES (1000,'a',1),(2000,'b',2),(3000,'a',3);
SELECT h AS hostname, sum(v) AS total FROM gap GROUP BY hostname ORDER BY hostname;
SELECT h, sum(v) FROM gap GROUP BY 1 ORDER BY 1;
SELECT h, sum(v) AS total FROM gap GROUP BY h ORDER BY total DESC;
