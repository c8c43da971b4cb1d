CREATE TABLE tb (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO tb (h, ts, v) VALUES ('a', 0, 1.0), ('a', 30000, 3.0), ('a', 60000, 5.0), ('b', 61000, 7.0);
SELECT date_trunc('minute', ts) AS m, max(v) FROM tb GROUP BY m ORDER BY m;
SELECT date_bin(INTERVAL '30 seconds', ts) AS b, count(*) FROM tb GROUP BY b ORDER BY b;
SELECT date_trunc('minute', ts) AS m, h, avg(v) FROM tb GROUP BY m, h ORDER BY m, h;
