CREATE TABLE tsq (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tsq VALUES (0,'a',0),(15000,'a',5),(30000,'a',20),(45000,'a',22),(60000,'a',50);
TQL EVAL (60, 60, '30s') max_over_time(sum(tsq)[1m:15s]);
TQL EVAL (60, 60, '30s') avg_over_time(rate(tsq[30s])[30s:15s]);
