CREATE TABLE tg (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tg VALUES (30000,'a',1),(30000,'b',0.5);
TQL EVAL (30, 30, '30s') sin(tg * 0) + cos(tg * 0);
TQL EVAL (30, 30, '30s') round(deg(tg), 0.001);
TQL EVAL (30, 30, '30s') round(atan(tg), 0.001);
TQL EVAL (30, 30, '30s') tg * 0 + pi();
TQL EVAL (30, 30, '30s') sort_by_label_desc(tg, 'h');
