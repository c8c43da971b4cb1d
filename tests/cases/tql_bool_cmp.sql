CREATE TABLE tbc (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tbc VALUES (30000,'a',1),(30000,'b',5),(30000,'c',9);
TQL EVAL (30, 30, '30s') tbc > 4;
TQL EVAL (30, 30, '30s') tbc > bool 4;
TQL EVAL (30, 30, '30s') tbc == 5;
TQL EVAL (30, 30, '30s') tbc != bool 5;
