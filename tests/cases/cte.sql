CREATE TABLE c1 (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO c1 (h, ts, v) VALUES ('a',1,1.0),('a',2,2.0),('b',1,3.0),('b',2,4.0);
WITH mx AS (SELECT h, max(v) AS mv FROM c1 GROUP BY h) SELECT h, mv FROM mx ORDER BY h;
WITH mx AS (SELECT h, max(v) AS mv FROM c1 GROUP BY h), big AS (SELECT h FROM mx WHERE mv > 2) SELECT count(*) AS c FROM big;
WITH x AS (SELECT 1 AS one) SELECT one + 1 AS two FROM x
