CREATE TABLE tm (h STRING, ts TIMESTAMP TIME INDEX, val DOUBLE, PRIMARY KEY (h));
INSERT INTO tm (h, ts, val) VALUES ('a',0,1.0),('a',15000,2.0),('a',30000,3.0),('b',0,10.0),('b',15000,20.0),('b',30000,30.0);
TQL EVAL (0, 30, '15s') tm;
TQL EVAL (30, 30, '15s') sum(tm);
TQL EVAL (30, 30, '15s') avg by (h) (tm)
