CREATE TABLE tsk (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tsk VALUES (30000,'a',5),(30000,'b',1),(30000,'c',9),(30000,'d',3);
TQL EVAL (30, 30, '30s') sort(tsk);
TQL EVAL (30, 30, '30s') sort_desc(tsk);
TQL EVAL (30, 30, '30s') topk(2, tsk);
TQL EVAL (30, 30, '30s') bottomk(2, tsk);
TQL EVAL (30, 30, '30s') quantile(0.5, tsk);
TQL EVAL (30, 30, '30s') count_values('band', tsk);
