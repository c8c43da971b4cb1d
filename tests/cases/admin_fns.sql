CREATE TABLE ad (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO ad (h, ts, v) VALUES ('a',1,1.0),('b',2,2.0);
ADMIN flush_table('ad');
SELECT count(*) FROM ad;
ADMIN compact_table('ad');
ADMIN gc(0);
SELECT count(*) FROM ad
