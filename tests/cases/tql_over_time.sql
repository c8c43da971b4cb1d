CREATE TABLE tot (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tot VALUES (0,'a',1),(15000,'a',3),(30000,'a',2),(45000,'a',8),(60000,'a',4);
TQL EVAL (60, 60, '30s') min_over_time(tot[1m]);
TQL EVAL (60, 60, '30s') max_over_time(tot[1m]);
TQL EVAL (60, 60, '30s') stddev_over_time(tot[1m]);
TQL EVAL (60, 60, '30s') quantile_over_time(0.5, tot[1m]);
TQL EVAL (60, 60, '30s') count_over_time(tot[1m]);
TQL EVAL (60, 60, '30s') last_over_time(tot[1m]);
