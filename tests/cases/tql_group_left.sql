CREATE TABLE tgl (ts TIMESTAMP TIME INDEX, job STRING, inst STRING, v DOUBLE, PRIMARY KEY (job, inst));
CREATE TABLE tglinfo (ts TIMESTAMP TIME INDEX, job STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tgl VALUES (30000,'api','i1',10),(30000,'api','i2',20),(30000,'db','i1',5);
INSERT INTO tglinfo VALUES (30000,'api',2),(30000,'db',4);
TQL EVAL (30, 30, '30s') tgl * on (job) group_left tglinfo;
TQL EVAL (30, 30, '30s') tgl / ignoring (inst) group_left tglinfo;
