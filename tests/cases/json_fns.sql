CREATE TABLE jf (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, doc STRING);
INSERT INTO jf VALUES (1000,'a','{"user":"u1","n":3,"ok":true}'),(2000,'b','{"user":"u2","n":7,"nested":{"x":1.5}}');
SELECT h, json_get_string(doc, '$.user') FROM jf ORDER BY h;
SELECT h, json_get_int(doc, '$.n') FROM jf ORDER BY h;
SELECT h, json_get_float(doc, '$.nested.x') FROM jf ORDER BY h;
SELECT h, json_path_exists(doc, '$.nested') FROM jf ORDER BY h;
