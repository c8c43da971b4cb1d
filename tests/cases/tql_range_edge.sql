CREATE TABLE tre (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tre VALUES (0,'a',10),(30000,'a',40);
TQL EVAL (0, 60, '30s') tre;
TQL EVAL (0, 60, '30s') rate(tre[30s]);
TQL EVAL (0, 60, '30s') increase(tre[1m]);
TQL EVAL (300, 330, '30s') tre;
