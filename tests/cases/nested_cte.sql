CREATE TABLE nc (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO nc VALUES (1000,'a',1),(2000,'b',2),(3000,'c',3),(4000,'a',4);
WITH s AS (SELECT h, sum(v) AS total FROM nc GROUP BY h), big AS (SELECT h, total FROM s WHERE total > 1) SELECT h, total FROM big ORDER BY total DESC;
WITH a AS (SELECT h, v FROM nc WHERE v > 1), b AS (SELECT h, v * 10 AS w FROM a) SELECT h, w FROM b ORDER BY w;
