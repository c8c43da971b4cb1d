CREATE TABLE he (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO he VALUES (1000,'a',1),(2000,'a',5),(3000,'b',2),(4000,'b',2),(5000,'c',9);
SELECT h, sum(v) FROM he GROUP BY h HAVING sum(v) > 4 ORDER BY h;
SELECT h, avg(v) FROM he GROUP BY h HAVING count(*) > 1 AND avg(v) < 4 ORDER BY h;
SELECT h, sum(v * 2) FROM he GROUP BY h HAVING sum(v * 2) >= 8 ORDER BY h;
