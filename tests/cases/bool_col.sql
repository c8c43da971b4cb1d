CREATE TABLE bc (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, ok BOOLEAN, v DOUBLE);
INSERT INTO bc VALUES (1000,'a',true,1),(2000,'b',false,2),(3000,'c',true,3);
SELECT h, ok FROM bc ORDER BY h;
SELECT count(*) FROM bc WHERE ok = true;
SELECT sum(v) FROM bc WHERE ok = false;
