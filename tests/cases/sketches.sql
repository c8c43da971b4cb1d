CREATE TABLE sk (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, u STRING, PRIMARY KEY (h)) WITH ('append_mode'='true');
INSERT INTO sk (h, ts, v, u) VALUES ('a',1,1.5,'u1'),('a',2,2.5,'u2'),('a',3,3.5,'u1'),('b',4,10.0,'u3'),('b',5,20.0,'u4');
SELECT approx_percentile(v, 0.5) AS med FROM sk;
SELECT h, approx_percentile(v, 1.0) AS mx FROM sk GROUP BY h ORDER BY h;
SELECT median(v) AS m FROM sk WHERE h = 'b';
SELECT hll_count(hll(u)) AS du FROM sk
