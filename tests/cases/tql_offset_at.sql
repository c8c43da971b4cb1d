CREATE TABLE toa (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO toa VALUES (0,'a',1),(15000,'a',2),(30000,'a',3),(45000,'a',4);
TQL EVAL (45, 45, '15s') toa;
TQL EVAL (45, 45, '15s') toa offset 15s;
TQL EVAL (45, 45, '15s') toa @ 15;
TQL EVAL (45, 45, '15s') toa offset 15s + toa;
