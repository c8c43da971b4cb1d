CREATE TABLE wf (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h));
INSERT INTO wf (h, ts, v) VALUES ('a',1,1.0),('a',2,2.0),('a',3,3.0),('b',1,10.0),('b',2,20.0);
SELECT h, ts, v, row_number() OVER (PARTITION BY h ORDER BY ts) AS rn FROM wf ORDER BY h, ts;
SELECT h, ts, sum(v) OVER (PARTITION BY h ORDER BY ts) AS run FROM wf ORDER BY h, ts;
SELECT h, ts, lag(v) OVER (PARTITION BY h ORDER BY ts) AS prev FROM wf ORDER BY h, ts
