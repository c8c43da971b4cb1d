CREATE TABLE ie (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO ie (h, ts, v) VALUES ('dup', 100, 1.0);
INSERT INTO ie (h, ts, v) VALUES ('dup', 100, 2.0);
SELECT h, ts, v FROM ie;
INSERT INTO ie (h, ts) VALUES ('noval', 200);
SELECT h, ts, v FROM ie ORDER BY ts;
INSERT INTO ie (ts, v, h) VALUES (300, 3.0, 'reorder');
SELECT h, v FROM ie WHERE ts = 300
