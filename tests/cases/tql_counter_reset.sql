CREATE TABLE tcr (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tcr VALUES (0,'a',100),(15000,'a',200),(30000,'a',10),(45000,'a',110),(60000,'a',210);
TQL EVAL (60, 60, '60s') rate(tcr[1m]);
TQL EVAL (60, 60, '60s') increase(tcr[1m]);
TQL EVAL (60, 60, '60s') resets(tcr[1m]);
