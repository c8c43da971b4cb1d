CREATE TABLE tqa (h STRING, ts TIMESTAMP TIME INDEX, val DOUBLE, PRIMARY KEY (h));
INSERT INTO tqa (h, ts, val) VALUES ('x',0,1.0),('x',15000,3.0),('y',0,2.0),('y',15000,6.0);
TQL EVAL (15, 15, '15s') sum(tqa);
TQL EVAL (15, 15, '15s') max(tqa);
TQL EVAL (15, 15, '15s') count(tqa);
TQL EVAL (15, 15, '15s') topk(1, tqa)
