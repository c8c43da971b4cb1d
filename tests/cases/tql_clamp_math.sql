CREATE TABLE tcm (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tcm VALUES (30000,'a',-2),(30000,'b',4),(30000,'c',9);
TQL EVAL (30, 30, '30s') clamp(tcm, 0, 5);
TQL EVAL (30, 30, '30s') clamp_min(tcm, 0);
TQL EVAL (30, 30, '30s') clamp_max(tcm, 5);
TQL EVAL (30, 30, '30s') abs(tcm);
TQL EVAL (30, 30, '30s') sgn(tcm);
TQL EVAL (30, 30, '30s') sqrt(clamp_min(tcm, 0));
TQL EVAL (30, 30, '30s') round(tcm / 4, 0.5);
