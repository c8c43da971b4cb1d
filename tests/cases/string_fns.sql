CREATE TABLE sf (h STRING, ts TIMESTAMP TIME INDEX, msg STRING, PRIMARY KEY (h));
INSERT INTO sf (h, ts, msg) VALUES ('a', 1, 'Hello World'), ('b', 2, 'greptime DB');
SELECT h, upper(msg) AS u FROM sf ORDER BY h;
SELECT h, lower(msg) AS l FROM sf ORDER BY h;
SELECT h, length(msg) AS n FROM sf ORDER BY h;
SELECT h, substr(msg, 1, 5) AS s FROM sf ORDER BY h;
SELECT h, concat(h, '-', msg) AS c FROM sf ORDER BY h;
SELECT h, replace(msg, 'o', '0') AS r FROM sf ORDER BY h;
SELECT h, trim('  pad  ') AS t FROM sf WHERE h = 'a'
