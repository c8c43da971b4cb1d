CREATE TABLE sm (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, s STRING);
INSERT INTO sm VALUES (1000,'a','Hello World'),(2000,'b','  pad  ');
SELECT h, upper(s), lower(s) FROM sm ORDER BY h;
SELECT h, length(s), char_length(s) FROM sm ORDER BY h;
SELECT h, substr(s, 1, 5) FROM sm ORDER BY h;
SELECT h, replace(s, 'l', 'L') FROM sm ORDER BY h;
SELECT h, concat(s, '!') FROM sm ORDER BY h;
SELECT h, trim(s) FROM sm ORDER BY h;
SELECT h, nullif(s, 'Hello World') FROM sm ORDER BY h;
