CREATE TABLE tab (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tab VALUES (0,'a',1),(30000,'a',2);
TQL EVAL (30, 30, '30s') absent(tab{h='zzz'});
TQL EVAL (30, 30, '30s') absent(tab{h='a'});
TQL EVAL (30, 30, '30s') absent_over_time(tab{h='zzz'}[1m]);
TQL EVAL (30, 30, '30s') absent_over_time(tab{h='a'}[1m]);
