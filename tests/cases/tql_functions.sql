CREATE TABLE tf (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tf VALUES (0,'a',0),(15000,'a',10),(30000,'a',20),(45000,'a',25),(60000,'a',50),(0,'b',5),(15000,'b',5),(30000,'b',2),(45000,'b',8),(60000,'b',8);
TQL EVAL (60, 60, '30s') deriv(tf[1m]);
TQL EVAL (60, 60, '30s') changes(tf[1m]);
TQL EVAL (60, 60, '30s') resets(tf[1m]);
TQL EVAL (60, 60, '30s') idelta(tf[1m]);
TQL EVAL (60, 60, '30s') predict_linear(tf[1m], 30);
