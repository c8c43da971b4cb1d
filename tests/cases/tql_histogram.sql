CREATE TABLE hq (ts TIMESTAMP TIME INDEX, le STRING, job STRING, v DOUBLE, PRIMARY KEY (job, le));
INSERT INTO hq VALUES (30000,'0.1','api',10),(30000,'0.5','api',30),(30000,'1','api',45),(30000,'+Inf','api',50);
INSERT INTO hq VALUES (30000,'0.1','db',5),(30000,'0.5','db',5),(30000,'1','db',20),(30000,'+Inf','db',20);
TQL EVAL (30, 30, '30s') histogram_quantile(0.5, hq);
TQL EVAL (30, 30, '30s') histogram_quantile(0.9, hq);
TQL EVAL (30, 30, '30s') histogram_quantile(0.99, sum by (le) (hq));
