CREATE TABLE rb (h STRING, r STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h, r));
INSERT INTO rb (h, r, ts, v) VALUES ('a','e',0,1.0),('a','e',10000,2.0),('b','w',0,5.0),('b','w',10000,6.0);
SELECT ts, h, sum(v) RANGE '10s' AS s FROM rb ALIGN '10s' BY (h) ORDER BY h, ts;
SELECT ts, max(v) RANGE '20s' AS m FROM rb ALIGN '10s' BY () ORDER BY ts
