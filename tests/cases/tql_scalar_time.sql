CREATE TABLE tst (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tst VALUES (0,'a',1),(15000,'a',3),(30000,'a',-2);
TQL EVAL (0, 30, '15s') time();
TQL EVAL (30, 30, '30s') tst * 0 + time();
TQL EVAL (30, 30, '30s') scalar(tst{h='a'}) * 2;
TQL EVAL (30, 30, '30s') vector(42);
TQL EVAL (30, 30, '30s') timestamp(tst) - time();
