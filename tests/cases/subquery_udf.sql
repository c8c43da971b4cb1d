CREATE TABLE su (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, doc JSON, PRIMARY KEY (h));
INSERT INTO su (h, ts, v, doc) VALUES ('a', 1000, 1.0, '{"k": 1}'), ('b', 2000, 5.0, '{"k": 2}'), ('c', 3000, 9.0, '{"nested": {"x": "y"}}');
SELECT h FROM su WHERE v > (SELECT avg(v) FROM su) ORDER BY h;
SELECT h, v FROM su WHERE h IN (SELECT h FROM su WHERE v >= 5.0) ORDER BY h;
SELECT h, v - (SELECT min(v) FROM su) AS d FROM su ORDER BY h;
SELECT upper(h) AS u, length(h) AS l FROM su ORDER BY h;
SELECT json_get_int(doc, 'k') AS k FROM su ORDER BY ts;
SELECT json_get_string(doc, 'nested.x') AS nx FROM su WHERE h = 'c';
SHOW DATABASES;
