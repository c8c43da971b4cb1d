CREATE TABLE dmr (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO dmr (h, ts, v) VALUES ('a',1,1.0),('a',2,2.0),('b',1,3.0);
DELETE FROM dmr WHERE h = 'a' AND ts = 1;
SELECT h, ts, v FROM dmr ORDER BY h, ts;
DELETE FROM dmr;
SELECT count(*) FROM dmr
