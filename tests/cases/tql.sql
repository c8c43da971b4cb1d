CREATE TABLE m1 (job STRING, ts TIMESTAMP TIME INDEX, greptime_value DOUBLE, PRIMARY KEY (job));
INSERT INTO m1 (job, ts, greptime_value) VALUES ('a', 0, 0.0), ('a', 30000, 30.0), ('a', 60000, 60.0), ('b', 60000, 5.0);
TQL EVAL (60, 60, '30s') m1;
TQL EVAL (60, 60, '30s') sum(m1);
TQL EVAL (60, 60, '30s') rate(m1{job="a"}[1m]);
